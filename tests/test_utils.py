"""Misc utils: memory probe string, Chinese char tokenization."""
from fengshen_amd.utils.utils import chinese_char_tokenize, report_memory


def test_report_memory_cpu_safe():
    s = report_memory("probe")
    assert "probe" in s and "memory" in s


def test_chinese_char_tokenize():
    out = chinese_char_tokenize("abc中文def")
    # CJK chars get space-separated; latin runs preserved
    assert "中" in out.split() and "文" in out.split()
    assert "abc" in out
    # idempotent-ish: already separated text keeps tokens
    again = chinese_char_tokenize(out)
    assert "中" in again.split()

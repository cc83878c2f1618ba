"""CLUE-1.1 task data -> UniMC unified format converters
(reference examples/clue1.1/data_preprocessing/*_preprocessing.py).

Each converter maps a CLUE jsonl line to
{"texta", "textb", "question", "choice", "answer", "label", "id"} —
the UniMC label-as-option input (clue1.1/solution/clue_unimc.py recipe).

Usage: python clue2unidata.py --task tnews --input train.json \
       --output tnews_train.jsonl
"""
from __future__ import annotations

import argparse
import json

TNEWS_LABELS = {
    "news_story": "故事", "news_culture": "文化",
    "news_entertainment": "娱乐", "news_sports": "体育",
    "news_finance": "财经", "news_house": "房产", "news_car": "汽车",
    "news_edu": "教育", "news_tech": "科技", "news_military": "军事",
    "news_travel": "旅游", "news_world": "国际", "news_stock": "股票",
    "news_agriculture": "农业", "news_game": "电竞",
}

IFLYTEK_QUESTION = "下面软件属于哪一个类别？"


def convert_tnews(data: dict) -> dict:
    choice = list(TNEWS_LABELS.values())
    answer = TNEWS_LABELS.get(data.get("label_desc", ""), "")
    return {
        "texta": data["sentence"], "textb": "",
        "question": "下面新闻属于哪一个类别？", "choice": choice,
        "answer": answer,
        "label": choice.index(answer) if answer else 0,
        "id": data.get("id", 0),
    }


def convert_afqmc(data: dict) -> dict:
    choice = ["不同", "相同"]
    label = int(data["label"]) if "label" in data else 0
    return {
        "texta": data["sentence1"], "textb": data["sentence2"],
        "question": "下面两个句子的语义是相同还是不同？", "choice": choice,
        "answer": choice[label] if "label" in data else "",
        "label": label, "id": data.get("id", 0),
    }


def convert_ocnli(data: dict) -> dict:
    mapping = {"entailment": "蕴含", "contradiction": "矛盾",
               "neutral": "中立"}
    choice = list(mapping.values())
    answer = mapping.get(data.get("label", ""), "")
    return {
        "texta": data["sentence1"], "textb": data["sentence2"],
        "question": "前提和假设是什么关系？", "choice": choice,
        "answer": answer,
        "label": choice.index(answer) if answer else 0,
        "id": data.get("id", 0),
    }


convert_cmnli = convert_ocnli


def convert_iflytek(data: dict) -> dict:
    # iflytek carries its label inventory in the data: use label_des
    choice = data.get("_choices") or []
    answer = data.get("label_des", "")
    return {
        "texta": data["sentence"], "textb": "",
        "question": IFLYTEK_QUESTION, "choice": choice,
        "answer": answer,
        "label": choice.index(answer) if answer in choice else 0,
        "id": data.get("id", 0),
    }


def convert_csl(data: dict) -> dict:
    choice = ["不是", "是"]
    label = int(data["label"]) if "label" in data else 0
    return {
        "texta": "；".join(data.get("keyword", [])),
        "textb": data["abst"],
        "question": "摘要的关键词是否全部为真实关键词？",
        "choice": choice,
        "answer": choice[label] if "label" in data else "",
        "label": label, "id": data.get("id", 0),
    }


def convert_wsc(data: dict) -> dict:
    target = data["target"]
    choice = ["指代", "不指代"]
    label = 0 if data.get("label", "true") == "true" else 1
    q = (f"句子中的代词“{target['span2_text']}”是否指代"
         f"“{target['span1_text']}”？")
    return {
        "texta": data["text"], "textb": "", "question": q,
        "choice": choice,
        "answer": choice[label] if "label" in data else "",
        "label": label, "id": data.get("id", 0),
    }


CONVERTERS = {
    "tnews": convert_tnews, "afqmc": convert_afqmc,
    "ocnli": convert_ocnli, "cmnli": convert_cmnli,
    "iflytek": convert_iflytek, "csl": convert_csl, "wsc": convert_wsc,
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--task", required=True, choices=sorted(CONVERTERS))
    p.add_argument("--input", required=True)
    p.add_argument("--output", required=True)
    args = p.parse_args()
    conv = CONVERTERS[args.task]
    with open(args.input, encoding="utf8") as f, \
            open(args.output, "w", encoding="utf8") as out:
        for line in f:
            if line.strip():
                out.write(json.dumps(conv(json.loads(line)),
                                     ensure_ascii=False) + "\n")


if __name__ == "__main__":
    main()

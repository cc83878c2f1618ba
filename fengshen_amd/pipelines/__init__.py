"""Task pipelines (reference fengshen/pipelines): each Pipeline wires a
model + tokenizer + collator + Trainer; `__call__` predicts, `.train()`
fine-tunes."""

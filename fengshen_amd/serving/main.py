"""FastAPI model server — one pipeline per process.

Behavioral parity: reference fengshen/API/main.py:12-74 (JSON config ->
pipeline import by type -> POST endpoint wrapping pipeline(input_text),
CORS, logging) + utils.py:14-128 (pydantic request/config).
Run: python -m fengshen_amd.serving.main config.json
"""
from __future__ import annotations

import json
import logging
import sys
from importlib import import_module
from typing import Any, Optional

from pydantic import BaseModel

logger = logging.getLogger("fengshen_amd.serving")


class RequestData(BaseModel):
    input_text: str
    extra: Optional[dict] = None


class GenerateRequest(BaseModel):
    input_text: str
    max_new_tokens: int = 64
    do_sample: bool = False
    top_k: int = 0
    top_p: float = 1.0
    temperature: float = 1.0


class APIConfig(BaseModel):
    pipeline_type: str = "text_classification"
    model: Optional[str] = None
    host: str = "127.0.0.1"
    port: int = 8000
    allow_origins: list = ["*"]
    pipeline_kwargs: dict = {}
    log_file: Optional[str] = None  # file+console logging (ref API/utils.py:131-155)

    @classmethod
    def from_json(cls, path: str) -> "APIConfig":
        with open(path) as f:
            return cls(**json.load(f))


def build_app(config: APIConfig, pipeline=None):
    from fastapi import FastAPI
    from fastapi.middleware.cors import CORSMiddleware

    if pipeline is None:
        mod = import_module(
            f"fengshen_amd.pipelines.{config.pipeline_type}")
        from transformers import AutoTokenizer
        tokenizer = AutoTokenizer.from_pretrained(config.model) \
            if config.model else None
        pipeline = mod.Pipeline(model=config.model, tokenizer=tokenizer,
                                **config.pipeline_kwargs)

    if config.log_file:
        fh = logging.FileHandler(config.log_file)
        fh.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)s %(message)s"))
        logger.addHandler(fh)
        logger.setLevel(logging.INFO)

    app = FastAPI(title=f"fengshen_amd {config.pipeline_type}")
    app.add_middleware(
        CORSMiddleware, allow_origins=config.allow_origins,
        allow_credentials=True, allow_methods=["*"], allow_headers=["*"])

    @app.get("/health")
    def health():
        return {"status": "ok", "pipeline": config.pipeline_type}

    @app.post("/predict")
    def predict(req: RequestData) -> Any:
        logger.info("request: %.80s", req.input_text)
        result = pipeline(req.input_text)
        return {"result": result}

    if hasattr(pipeline, "generate"):
        # generation endpoint (e.g. a model wrapped with the hipGraph
        # GraphedDecoder for launch-free decode; ref alt serving path
        # examples/mt5_summary/fastapi_mt5_summary.py:28-40)
        @app.post("/generate")
        def generate(req: GenerateRequest) -> Any:
            logger.info("generate: %.80s", req.input_text)
            kw = {}
            if req.do_sample:
                kw = dict(do_sample=True, top_k=req.top_k,
                          top_p=req.top_p, temperature=req.temperature)
            out = pipeline.generate(req.input_text,
                                    max_new_tokens=req.max_new_tokens, **kw)
            return {"result": out}

    return app


def main():
    cfg_path = sys.argv[1] if len(sys.argv) > 1 else None
    config = APIConfig.from_json(cfg_path) if cfg_path else APIConfig()
    app = build_app(config)
    import uvicorn
    uvicorn.run(app, host=config.host, port=config.port)


if __name__ == "__main__":
    main()

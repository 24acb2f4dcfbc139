"""RAG eval-suite driver.

Reference parity: distllm/rag/evaluate.py:18-99
(RetrievalAugmentedGenerationConfig, EvalSuiteConfig, run_eval_suite).
"""

from __future__ import annotations

import argparse
import json
from pathlib import Path
from typing import Any

from pydantic import Field

from ..generate import get_generator
from ..utils import BaseConfig
from .response_synthesizer import RagGenerator
from .search import RetrieverConfig
from .tasks import get_task


class RetrievalAugmentedGenerationConfig(BaseConfig):
    generator_config: dict[str, Any]
    retriever_config: RetrieverConfig | None = None

    def get_rag_model(self) -> RagGenerator:
        generator = get_generator(dict(self.generator_config))
        retriever = (
            self.retriever_config.get_retriever()
            if self.retriever_config is not None
            else None
        )
        return RagGenerator(generator=generator, retriever=retriever)


class EvalSuiteConfig(BaseConfig):
    rag_configs: list[RetrievalAugmentedGenerationConfig]
    tasks: list[str] = Field(description="task names (see rag.tasks.TASKS)")
    download_dir: Path = Path("eval_data")
    output_path: Path = Path("eval_results.json")


def run_eval_suite(config: EvalSuiteConfig) -> list[dict[str, Any]]:
    results = []
    for rag_config in config.rag_configs:
        model = rag_config.get_rag_model()
        for task_name in config.tasks:
            task = get_task(task_name, config.download_dir)
            scores = task.evaluate(model)
            row = {
                "task": task_name,
                "generator": rag_config.generator_config.get("name"),
                "rag": rag_config.retriever_config is not None,
                **scores,
            }
            print(row)
            results.append(row)
    config.output_path.parent.mkdir(parents=True, exist_ok=True)
    config.output_path.write_text(json.dumps(results, indent=2))
    return results


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--config", type=Path, required=True)
    args = parser.parse_args()
    run_eval_suite(EvalSuiteConfig.from_yaml(args.config))

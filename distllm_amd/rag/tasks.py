"""RAG evaluation tasks (QA benchmarks).

Reference parity: distllm/rag/tasks/* (litqa, sciq, pubmedqa,
protein_function_qa, protein_interaction_qa). Same data formats,
download URLs and MC-question construction; download uses curl (this
environment is offline — pre-place the files under download_dir to run).

Deviation noted: the reference's compute_precision zips ground truths
against the *filtered* predictions (misaligned pairs,
rag/tasks/base.py:107-130); here abstained pairs are dropped together.
"""

from __future__ import annotations

import json
import random
import subprocess
from abc import ABC, abstractmethod
from pathlib import Path

from ..generate.prompts import get_prompt_template

ABSTAIN = "I cannot answer."


def curl_download(out_path: Path, url: str) -> None:
    if Path(out_path).exists():
        return
    subprocess.run(["curl", "-L", "-o", str(out_path), url], check=True)


class QuestionAnswerTask(ABC):
    task_name = ""
    download_url = ""

    def __init__(self, download_dir: Path):
        if not self.task_name:
            raise NotImplementedError("task_name must be set")
        self.prompt_template = get_prompt_template(
            {"name": "question_answer"}
        )
        self.download_dir = Path(download_dir) / self.task_name
        self.download_dir.mkdir(parents=True, exist_ok=True)
        self.data_file = self.download_dir / Path(self.download_url).name

    def download(self) -> None:
        curl_download(self.data_file, self.download_url)

    @abstractmethod
    def load_data(self) -> tuple[list[str], list[str]]:
        ...

    @staticmethod
    def _mc(question: str, options: list[str]) -> str:
        mark = "" if question.endswith("?") else "?"
        lines = [f"{question}{mark}", "Options:"]
        lines += [f"{i + 1}. {o}" for i, o in enumerate(options)]
        return "\n".join(lines) + "\n"

    def compute_accuracy(self, ground_truths, preds) -> float:
        if not ground_truths:
            return 0.0
        correct = sum(g == a for g, a in zip(ground_truths, preds))
        return correct / len(ground_truths)

    def compute_precision(self, ground_truths, preds) -> float:
        pairs = [(g, a) for g, a in zip(ground_truths, preds)
                 if a != ABSTAIN]
        if not pairs:
            return 0.0
        return sum(g == a for g, a in pairs) / len(pairs)

    def evaluate(self, generator) -> dict[str, float]:
        self.download()
        questions, ground_truths = self.load_data()
        preds = generator.generate(questions, self.prompt_template)
        return {
            "accuracy": self.compute_accuracy(ground_truths, preds),
            "precision": self.compute_precision(ground_truths, preds),
        }


class LitQATask(QuestionAnswerTask):
    task_name = "litqa"
    download_url = ("https://raw.githubusercontent.com/Future-House/"
                    "LitQA/main/litqa-v0.jsonl")

    def load_data(self):
        questions, answers = [], []
        for line in self.data_file.read_text().splitlines():
            if not line.strip():
                continue
            e = json.loads(line)
            ideal = e["ideal"].lower()
            distractors = [d.lower() for d in e["distractors"]]
            chosen = random.sample(distractors, min(3, len(distractors)))
            chosen += [""] * (3 - len(chosen))
            options = [ideal, *chosen]
            random.shuffle(options)
            questions.append(self._mc(e["question"], options))
            answers.append(ideal)
        return questions, answers


class SciQTask(QuestionAnswerTask):
    task_name = "sciq"
    download_url = ("https://raw.githubusercontent.com/ogkdmr/"
                    "sciqa_questions/main/test.json")

    def load_data(self):
        data = json.loads(self.data_file.read_text())
        questions, answers = [], []
        for e in data:
            options = [e["correct_answer"], e["distractor1"],
                       e["distractor2"], e["distractor3"]]
            questions.append(self._mc(e["question"], options))
            answers.append(e["correct_answer"].lower())
        return questions, answers


class PubmedQATask(QuestionAnswerTask):
    task_name = "pubmedqa"
    download_url = ("https://raw.githubusercontent.com/pubmedqa/"
                    "pubmedqa/master/data/ori_pqal.json")

    def load_data(self):
        data = json.loads(self.data_file.read_text())
        questions, answers = [], []
        for _k, e in data.items():
            q = e["QUESTION"]
            mark = "" if q.endswith("?") else "?"
            ctx = "\n".join(e["CONTEXTS"])
            mc = (f"Most relevant context:\n{ctx}\n{q}{mark}\n"
                  "Options:\n1. yes\n2. no\n3. maybe\n")
            questions.append(mc)
            answers.append(e["final_decision"].lower())
        return questions, answers


class ProteinFunctionQATask(QuestionAnswerTask):
    task_name = "protein_function_qa"
    download_url = ("https://raw.githubusercontent.com/ramanathanlab/"
                    "AmpQA/main/FunctionQA.jsonl")
    max_question_words = 200

    def load_data(self):
        questions, answers = [], []
        for line in self.data_file.read_text().splitlines():
            if not line.strip():
                continue
            e = json.loads(line)
            q = e.get("question", "")
            if len(q.split()) > self.max_question_words:
                continue
            options = [str(o) for o in e.get("options", [])]
            if options:
                questions.append(self._mc(q, options))
            else:
                questions.append(q)
            answers.append(str(e.get("answer", "")).lower())
        return questions, answers


class ProteinInteractionQATask(QuestionAnswerTask):
    task_name = "protein_interaction_qa"
    download_url = ("https://raw.githubusercontent.com/ramanathanlab/"
                    "AmpQA/main/interactionQA.json")

    def load_data(self):
        data = json.loads(self.data_file.read_text())
        entries = data if isinstance(data, list) else [
            e for v in data.values() for e in (v if isinstance(v, list)
                                               else [v])
        ]
        questions, answers = [], []
        for e in entries:
            q = e.get("question", "")
            options = [str(o) for o in e.get("options", [])]
            questions.append(self._mc(q, options) if options else q)
            answers.append(str(e.get("answer", "")).lower())
        return questions, answers


TASKS: dict[str, type[QuestionAnswerTask]] = {
    "litqa": LitQATask,
    "sciq": SciQTask,
    "pubmedqa": PubmedQATask,
    "protein_function_qa": ProteinFunctionQATask,
    "protein_interaction_qa": ProteinInteractionQATask,
}


def get_task(name: str, download_dir: Path) -> QuestionAnswerTask:
    if name not in TASKS:
        raise ValueError(f"unknown task {name!r}; options: {sorted(TASKS)}")
    return TASKS[name](download_dir)

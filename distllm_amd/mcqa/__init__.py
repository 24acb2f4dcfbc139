from .harness import MCQAConfig, run_mcqa  # noqa: F401

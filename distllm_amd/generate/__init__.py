"""Generate subsystem: readers, prompts, generators, writers."""

from .generators import GeneratorConfigs, get_generator  # noqa: F401
from .prompts import PromptConfigs, get_prompt_template  # noqa: F401
from .readers import ReaderConfigs, get_reader  # noqa: F401
from .writers import GenWriterConfigs, get_generate_writer  # noqa: F401

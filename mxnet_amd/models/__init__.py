"""Model families beyond the vision model zoo (BERT, language models)."""
from . import bert  # noqa: F401

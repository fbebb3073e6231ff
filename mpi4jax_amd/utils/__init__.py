from .tokens import NOTSET, Token, raise_if_token_is_set  # noqa: F401
from .validation import enforce_types  # noqa: F401
from .status import Status, ANY_SOURCE, ANY_TAG  # noqa: F401

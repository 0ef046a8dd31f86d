from .token import CachedToken, TokenError  # noqa: F401
from .cm import FTICMClient  # noqa: F401
from .fm import FTIFMClient  # noqa: F401

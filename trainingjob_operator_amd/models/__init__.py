
# importing the package registers every model family in config.CONFIGS
from . import moe_llama  # noqa: E402,F401

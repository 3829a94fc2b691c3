from .containers import AttrDict
from .seed import fix_seed
from . import rotate

__all__ = ["AttrDict", "fix_seed", "rotate"]

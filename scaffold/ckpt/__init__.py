from .layout import latest_checkpoint, read_manifest, LAYOUT_VERSION
from .sharded import save_sharded, load_sharded

__all__ = ["latest_checkpoint", "read_manifest", "LAYOUT_VERSION", "save_sharded", "load_sharded"]

from .meta_allocator import (init_meta_allocator, load_allocator_ext,  # noqa: F401
                             allocator_installed)

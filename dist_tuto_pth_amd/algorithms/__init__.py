from .ring import ring_all_reduce, chunked_ring_all_reduce, gather_to_root  # noqa: F401

from .collectives import gather_embeddings, gather_ids, sync_sparse_grads, world_info

__all__ = ["gather_embeddings", "gather_ids", "sync_sparse_grads", "world_info"]

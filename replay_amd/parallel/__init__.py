from .collectives import gather_embeddings, gather_ids, world_info

__all__ = ["gather_embeddings", "gather_ids", "world_info"]

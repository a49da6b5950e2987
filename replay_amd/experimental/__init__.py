"""Experimental namespace (reference replay/experimental — layer 12)."""

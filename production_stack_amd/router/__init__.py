"""Router package (filled in router milestone)."""

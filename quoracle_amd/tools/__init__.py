"""Dev tooling (the reference's mix tasks)."""

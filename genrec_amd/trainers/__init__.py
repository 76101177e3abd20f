"""Trainer entry points.

Each trainer module registers its own gin-configurable ``train`` — import
only the trainer you run (as the reference does via per-script configs);
importing several in one process re-registers the name, last one wins.
"""

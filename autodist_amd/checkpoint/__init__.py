"""Checkpointing (reference: autodist/checkpoint/)."""
from autodist_amd.checkpoint.saved_model_builder import SavedModelBuilder
from autodist_amd.checkpoint.saver import Saver

__all__ = ["Saver", "SavedModelBuilder"]

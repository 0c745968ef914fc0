"""Workflow runtime: train / deploy / batch-predict / eval executables."""

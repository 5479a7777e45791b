"""Default export generator (reference default_export_generator.py:33-133)."""

from __future__ import annotations

from tensor2robot_amd import gin
from tensor2robot_amd.export_generators import abstract_export_generator


@gin.configurable
class DefaultExportGenerator(abstract_export_generator.AbstractExportGenerator):
  """Numpy + tf_example receiver export (the standard servable)."""

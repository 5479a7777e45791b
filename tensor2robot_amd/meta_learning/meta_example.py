"""MetaExample packing: merge per-episode examples into one record.

Reference `meta_learning/meta_example.py:27-65`: make_meta_example packs
condition/inference episode tf.Examples into a single Example (or
SequenceExample) whose keys carry 'condition_ep{i}/' / 'inference_ep{i}/'
prefixes; the FixedLenMetaExamplePreprocessor's exploded spec
(preprocessors.create_metaexample_spec) parses them back.

Examples here are the native dict representation used by
tensor2robot_amd.data.example (feature dict, or (context, feature_lists)
tuple for sequence examples).
"""

from __future__ import annotations

from typing import Dict, List, Sequence, Tuple, Union

ExampleDict = Dict[str, object]
SequenceExampleTuple = Tuple[ExampleDict, Dict[str, Sequence[object]]]


def append_example(meta: ExampleDict, ep_example: ExampleDict,
                   prefix: str) -> None:
  """Adds an episode Example's features under prefix/ (reference :46-52)."""
  for key, value in ep_example.items():
    meta[f"{prefix}/{key}"] = value


def append_sequence_example(meta: SequenceExampleTuple,
                            ep_example: SequenceExampleTuple,
                            prefix: str) -> None:
  """Adds a SequenceExample's context + feature lists (reference :54-65)."""
  context, feature_lists = meta
  ep_context, ep_lists = ep_example
  for key, value in ep_context.items():
    context[f"{prefix}/{key}"] = value
  for key, value in ep_lists.items():
    feature_lists[f"{prefix}/{key}"] = value


def make_meta_example(
    condition_examples: List[Union[ExampleDict, SequenceExampleTuple]],
    inference_examples: List[Union[ExampleDict, SequenceExampleTuple]]):
  """Creates a single MetaExample (reference :27-43)."""
  first = condition_examples[0]
  is_sequence = isinstance(first, tuple)
  if is_sequence:
    meta: SequenceExampleTuple = ({}, {})
    append_fn = append_sequence_example
  else:
    meta = {}
    append_fn = append_example
  for i, ex in enumerate(condition_examples):
    append_fn(meta, ex, f"condition_ep{i}")
  for i, ex in enumerate(inference_examples):
    append_fn(meta, ex, f"inference_ep{i}")
  return meta

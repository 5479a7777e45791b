"""Run-mode constants (the Estimator ModeKeys analog)."""

TRAIN = "train"
EVAL = "eval"
PREDICT = "predict"

ALL_MODES = (TRAIN, EVAL, PREDICT)


def validate(mode):
  if mode not in ALL_MODES:
    raise ValueError(f"Unknown mode {mode!r}; expected one of {ALL_MODES}")
  return mode

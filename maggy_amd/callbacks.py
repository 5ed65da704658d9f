"""Training-loop callbacks that report metrics to the driver.

Parity: /root/reference/maggy/callbacks.py:20-66 — the reference ships
Keras ``on_batch_end``/``on_epoch_end`` callbacks; this framework is
PyTorch-only, so the same capability is a pair of small helpers for plain
torch training loops, plus duck-typed Keras-style classes for users porting
reference code.
"""


class BatchEnd:
    """Report a chosen metric every batch: ``cb = BatchEnd(reporter,
    "loss"); cb(step, {"loss": ...})``."""

    def __init__(self, reporter, metric="loss"):
        self.reporter = reporter
        self.metric = metric

    def __call__(self, step, logs):
        if logs and self.metric in logs:
            self.reporter.broadcast(logs[self.metric], step)


class EpochEnd:
    """Report a chosen metric once per epoch."""

    def __init__(self, reporter, metric="val_loss"):
        self.reporter = reporter
        self.metric = metric

    def __call__(self, epoch, logs):
        if logs and self.metric in logs:
            self.reporter.broadcast(logs[self.metric], epoch)


# Keras-style duck-typed names kept for reference-code ports
class KerasBatchEnd(BatchEnd):
    def on_batch_end(self, batch, logs=None):
        self(batch, logs or {})


class KerasEpochEnd(EpochEnd):
    def on_epoch_end(self, epoch, logs=None):
        self(epoch, logs or {})

"""Trainer callbacks.

Reference parity: pipegoose/trainer/callback.py:9-13 declared two events and
no dispatch; here the full event set is real and the Trainer fires them.
Callbacks are ordered by ``order`` (same convention as the pipeline job
callbacks, reference _job/callback.py:5-30).
"""


class Callback:
    order: int = 0

    def on_fit_start(self, trainer):        ...
    def on_fit_end(self, trainer):          ...
    def on_epoch_start(self, trainer):      ...
    def on_epoch_end(self, trainer):        ...
    def on_step_start(self, trainer):       ...
    def on_step_end(self, trainer, loss):   ...
    def on_eval_start(self, trainer):       ...
    def on_eval_end(self, trainer, metrics): ...


class CallbackList:
    def __init__(self, callbacks):
        self.callbacks = sorted(callbacks or [], key=lambda c: c.order)

    def fire(self, event: str, *args):
        for cb in self.callbacks:
            getattr(cb, event)(*args)

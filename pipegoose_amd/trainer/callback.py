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


class CheckpointCallback(Callback):
    """Periodic checkpoint: weights (async D2H+write) + optimizer/scheduler/
    step state in the reference's shard layout every ``every_steps`` steps
    and at fit end (nn/utils.py)."""

    def __init__(self, ckpt_dir: str, every_steps: int = 0):
        self.ckpt_dir = ckpt_dir
        self.every_steps = every_steps

    def _save(self, trainer):
        from pipegoose_amd.nn.utils import save_pretrained, save_training_state
        save_pretrained(trainer.model, self.ckpt_dir,
                        parallel_context=trainer.ctx, async_save=True)
        save_training_state(trainer.optimizer, self.ckpt_dir,
                            parallel_context=trainer.ctx,
                            step=trainer.state.global_step,
                            lr_scheduler=trainer.lr_scheduler)

    def on_step_end(self, trainer, loss):
        if self.every_steps and trainer.state.global_step % self.every_steps == 0:
            self._save(trainer)

    def on_fit_end(self, trainer):
        from pipegoose_amd.nn.utils import wait_for_async_saves
        self._save(trainer)
        wait_for_async_saves()

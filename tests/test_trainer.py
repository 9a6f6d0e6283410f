"""Trainer loop + callbacks + logger (real implementations of the reference's
trainer stubs, pipegoose/trainer/trainer.py:29-35)."""
import torch

from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
from pipegoose_amd.testing.utils import init_parallel_context, spawn
from pipegoose_amd.trainer import (Callback, DistributedLogger, ThroughputMeter,
                                   Trainer, TrainerStage)


class _Recorder(Callback):
    def __init__(self):
        self.events = []

    def on_fit_start(self, trainer):
        self.events.append("fit_start")

    def on_epoch_start(self, trainer):
        self.events.append("epoch_start")

    def on_step_end(self, trainer, loss):
        self.events.append(("step", trainer.state.global_step, loss))

    def on_fit_end(self, trainer):
        self.events.append("fit_end")


def _batches(n, batch=2, seq=16, vocab=256):
    torch.manual_seed(0)
    for _ in range(n):
        ids = torch.randint(0, vocab, (batch, seq))
        yield {"input_ids": ids, "labels": ids}


def _run_trainer_loop(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(5)
    model = BloomForCausalLM(bloom_tiny(), ctx)
    optim = torch.optim.SGD(model.parameters(), lr=0.05)
    rec = _Recorder()
    trainer = Trainer(model, optim, parallel_context=ctx, callbacks=[rec],
                      max_grad_norm=1.0, log_interval=0)
    state = trainer.fit(list(_batches(8)), epochs=1)

    assert state.stage == TrainerStage.FINISHED
    assert state.global_step == 8
    assert state.tokens_seen == 8 * 2 * 16
    assert rec.events[0] == "fit_start" and rec.events[-1] == "fit_end"
    steps = [e for e in rec.events if isinstance(e, tuple)]
    assert len(steps) == 8
    # training on repeated tiny data must reduce the loss
    assert steps[-1][2] < steps[0][2]

    metrics = trainer.evaluate(list(_batches(2)))
    assert "eval_loss" in metrics and metrics["eval_loss"] > 0
    ctx.destroy()


def test_trainer_fit_and_eval():
    spawn(_run_trainer_loop, world_size=1)


def test_throughput_meter():
    m = ThroughputMeter(window=4)
    for _ in range(6):
        m.update(100)
    assert m.tokens_per_sec > 0
    assert m.ms_per_step >= 0


def test_logger_rank_prefix(capsys):
    log = DistributedLogger("t_logger", parallel_context=None)
    log.info("hello")  # no context → always logs, no crash


def _run_trainer_with_pp2(rank, world_size, port):
    """Trainer driving a pipeline-parallel model (the examples path)."""
    from pipegoose_amd.models.bloom import (BloomForCausalLM, bloom_tiny,
                                            make_causal_lm_loss)
    from pipegoose_amd.nn.pipeline_parallel import PipelineParallel
    ctx = init_parallel_context(rank, world_size, port,
                                pipeline_parallel_size=2)
    torch.manual_seed(6)
    model = BloomForCausalLM(bloom_tiny(), ctx)
    model = PipelineParallel(model, ctx, n_microbatches=2,
                             loss_fn=make_causal_lm_loss(ctx)).parallelize()
    optim = torch.optim.SGD(model.parameters(), lr=0.01)
    trainer = Trainer(model, optim, parallel_context=ctx, log_interval=0)

    torch.manual_seed(7)
    batches = [{"input_ids": torch.randint(0, 256, (4, 16)),
                "labels": torch.randint(0, 256, (4, 16))} for _ in range(3)]
    # PP: loss returned only on the last stage; trainer must tolerate None
    state = trainer.fit(batches, epochs=1)
    assert state.global_step == 3
    ctx.destroy()


def test_trainer_with_pipeline_parallel():
    spawn(_run_trainer_with_pp2, world_size=2)

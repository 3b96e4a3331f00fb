"""Trainer metric publishing: the "<tag> last" loss must be the true
last-batch loss, not the window average again (reference trainer.py
publishes the last batch loss)."""

import torch

from modalities_amd.batch import DatasetBatch
from modalities_amd.logging_broker.broker import (MessageBroker, MessagePublisher,
                                                  MessageTypes)
from modalities_amd.loss_functions import CLMCrossEntropyLoss
from modalities_amd.training.trainer import Trainer


class _CaptureSubscriber:
    def __init__(self):
        self.messages = []

    def consume_message(self, message):
        self.messages.append(message)

    def consume_dict(self, message_dict):
        pass


class _TinyModel(torch.nn.Module):
    def __init__(self, vocab=17):
        super().__init__()
        self.emb = torch.nn.Embedding(vocab, 8)
        self.head = torch.nn.Linear(8, vocab)

    def forward(self, inputs):
        return {"logits": self.head(self.emb(inputs["input_ids"]))}


def _batches(n, vocab=17, seed=0):
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n):
        x = torch.randint(0, vocab, (2, 6), generator=g)
        out.append(DatasetBatch(samples={"input_ids": x[:, :-1]},
                                targets={"target_ids": x[:, 1:]}))
    return out


def test_last_loss_is_last_batch_not_window_average():
    torch.manual_seed(7)
    model = _TinyModel()
    opt = torch.optim.SGD(model.parameters(), lr=0.5)  # big lr -> losses move
    broker = MessageBroker()
    cap = _CaptureSubscriber()
    broker.add_subscriber(MessageTypes.EVALUATION_RESULT, cap)
    pub = MessagePublisher(broker, global_rank=0, local_rank=0)

    n_steps = 4
    trainer = Trainer(
        global_rank=0, progress_publisher=pub, evaluation_result_publisher=pub,
        gradient_acc_steps=1, global_num_tokens_per_train_step=10,
        num_seen_train_steps=0, global_num_seen_tokens=0,
        num_target_steps=n_steps, num_target_tokens=10 * n_steps,
        training_log_interval_in_steps=n_steps)  # one window over all steps

    loss_fun = CLMCrossEntropyLoss(target_key="target_ids",
                                   prediction_key="logits")
    trainer.train(model, _batches(n_steps), opt, None, loss_fun)

    evals = [m.payload for m in cap.messages
             if getattr(m.payload, "dataloader_tag", None) == "train"]
    assert evals, "no train metrics published"
    losses = evals[-1].losses
    avg = losses[f"{loss_fun.tag} average"].value
    last = losses[f"{loss_fun.tag} last"].value
    # with a large LR over 4 batches the last loss differs from the average
    assert abs(float(avg) - float(last)) > 1e-4, (avg, last)

    # and "last" matches an independent recompute of the final batch's loss
    torch.manual_seed(7)
    model2 = _TinyModel()
    opt2 = torch.optim.SGD(model2.parameters(), lr=0.5)
    seen = []
    for batch in _batches(n_steps):
        out = model2(batch.samples)
        loss = torch.nn.functional.cross_entropy(
            out["logits"].reshape(-1, 17), batch.targets["target_ids"].reshape(-1))
        seen.append(loss.item())
        loss.backward()
        opt2.step()
        opt2.zero_grad()
    assert float(last) == torch.tensor(seen[-1]).item()

"""Sequential (reference-faithful) local training engine: one client at a
time (reference: src/train_classifier_fed.py:106-121, 184-210).  This is the
semantics oracle; the MI355X fast path lives in fed/batched.py.
"""
import torch

from ..models import make_model
from ..utils import make_optimizer, collate, to_device
from ..metrics import Metric


class SequentialClientTrainer:
    """Trains one client's slice for cfg['num_epochs']['local'] epochs of
    momentum-SGD with global-norm grad clipping at 1.0."""

    def __init__(self, cfg):
        self.cfg = cfg
        self._model_cache = {}

    def _model_for(self, rate, device):
        key = (rate, str(device))
        if key not in self._model_cache:
            self._model_cache[key] = make_model(self.cfg, model_rate=rate).to(device)
        return self._model_cache[key]

    def train_clients(self, client_slots, user_idx, local_parameters,
                      model_rate, make_loader, label_split, lr, logger=None):
        """Train the given slots (indices into user_idx) one at a time.
        Returns [(slot, trained_state_dict), ...]."""
        out = []
        for m in client_slots:
            user = user_idx[m]
            loader = make_loader(user)
            trained = self.train_client(
                local_parameters[m], model_rate[user], loader,
                label_split[user], lr, logger)
            out.append((m, trained))
        return out

    def train_client(self, local_parameters, rate, data_loader, label_split,
                     lr, logger=None, metric=None, is_lm=False):
        cfg = self.cfg
        device = cfg['device']
        model = self._model_for(rate, device)
        model.load_state_dict(local_parameters)
        model.train(True)
        optimizer = make_optimizer(model, lr, cfg)
        metric = metric or Metric()
        for _ in range(cfg['num_epochs']['local']):
            for input in data_loader:
                input = collate(input)
                input_size = input['label'].size(0)
                input['label_split'] = torch.tensor(label_split)
                input = to_device(input, device)
                optimizer.zero_grad()
                output = model(input)
                output['loss'].backward()
                torch.nn.utils.clip_grad_norm_(model.parameters(), 1)
                optimizer.step()
                if logger is not None:
                    evaluation = metric.evaluate(
                        cfg['metric_name']['train']['Local'], input, output)
                    logger.append(evaluation, 'train', n=input_size)
        return {k: v.detach().clone() for k, v in model.state_dict().items()}

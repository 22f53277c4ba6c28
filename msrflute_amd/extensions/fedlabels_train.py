"""FedLabels semi-supervised client training.

Reference: core/trainer.py:503-619 (``run_train_epoch_sup``) plus the VAT
label-estimation helper (utils/utils.py:618-678).  Supervised warm-up on
the labeled shard, then (after the burn-out round) pseudo-label estimation
by comparing local vs. server logits and an unsupervised update of a
separate network trained with CE + KL-consistency + L2-to-initial losses.
"""

from __future__ import annotations

import copy
import random

import torch
from torch.utils.data import DataLoader, Subset

from ..utils import to_device


def get_label_VAT(local_logits, server_logits, thre, comp):
    """Estimate pseudo-labels by confidence/variance comparison
    (reference: utils/utils.py:618-678)."""
    bs = local_logits.shape[0]
    labels, idx, var = [], [], []
    server = local = 0
    ratio = 0

    if comp == "var":
        local_var = torch.var(local_logits, dim=1)
        server_var = torch.var(server_logits, dim=1)
        for i in range(bs):
            if local_var[i] >= server_var[i] and torch.max(local_logits[i]) > thre:
                labels.append(torch.argmax(local_logits[i]))
                idx.append(i)
                var.append(server_var[i] / local_var[i])
                local += 1
            if local_var[i] < server_var[i] and torch.max(server_logits[i]) > thre:
                labels.append(torch.argmax(server_logits[i]))
                idx.append(i)
                var.append(local_var[i] / server_var[i])
                server += 1
    elif comp == "ent":
        import scipy.stats as scipyst
        local_var = scipyst.entropy(local_logits.cpu(), axis=1) + 1e-5
        server_var = scipyst.entropy(server_logits.cpu(), axis=1) + 1e-5
        for i in range(bs):
            if 1 / local_var[i] >= 1 / server_var[i] and torch.max(local_logits[i]) > thre:
                labels.append(torch.argmax(local_logits[i]))
                idx.append(i)
                var.append(torch.tensor((1 / server_var[i]) / (1 / local_var[i])))
                local += 1
            if 1 / local_var[i] < 1 / server_var[i] and torch.max(server_logits[i]) > thre:
                labels.append(torch.argmax(server_logits[i]))
                idx.append(i)
                var.append(torch.tensor((1 / local_var[i]) / (1 / server_var[i])))
                server += 1

    if labels:
        labels = torch.stack(labels)
        var = torch.stack([torch.as_tensor(v) for v in var])
        ratio = server / (server + local)
    return labels, idx, var, ratio


def run_train_epoch_sup(trainer, desired_max_samples=None,
                        apply_privacy_metrics=False, algo_payload=None):
    """Returns (total_est_labels, train_loss, unsup_state_dict)."""
    sum_train_loss = 0.0
    num_samples = 0
    round_ = algo_payload["iter"]
    cfg = algo_payload["config"]
    trainer.reset_gradient_power()
    trainer.zero_grad()

    KL_pointLoss = torch.nn.KLDivLoss(reduction="none", log_target=True)
    MSELoss = torch.nn.MSELoss()
    LogSoftmax = torch.nn.LogSoftmax(dim=1)
    Softmax = torch.nn.Softmax(dim=1)
    loss_func = torch.nn.CrossEntropyLoss()
    initial_net = copy.deepcopy(trainer.model)

    normal_dataset, unsupdataset, unsupdataset_rand = algo_payload["data"]
    trainer.optimizer = torch.optim.SGD(trainer.model.parameters(), lr=0.003,
                                        momentum=0)

    # supervised warm-up (reference: trainer.py:535-546)
    loss = None
    for _ in range(int(cfg["train_ep"])):
        sup_train = DataLoader(normal_dataset, batch_size=64, shuffle=True)
        images, labels = next(iter(sup_train))
        trainer.zero_grad()
        labels = to_device(labels)
        log_probs = trainer.model(to_device(images))
        loss = loss_func(log_probs, labels)
        num_samples += len(labels)
        sum_train_loss += loss.item()
        loss.backward()
        trainer.optimizer.step()

    trainer.estimate_sufficient_stats()
    trainer.step += 1

    # unsupervised phase (reference: trainer.py:552-617)
    net = copy.deepcopy(initial_net)
    optimizer = torch.optim.SGD(net.parameters(), lr=cfg["eta"], momentum=0)
    total_est_labels = 0

    if round_ >= cfg["burnout_round"]:
        for _ in range(int(cfg["unsuptrain_ep"])):
            unl_bs = min(int(cfg["unl_bs"]), len(unsupdataset))
            if unl_bs == 0:
                break
            data_idx = random.sample(range(len(unsupdataset)), unl_bs)
            ldr = DataLoader(Subset(unsupdataset, indices=data_idx),
                             batch_size=cfg["bs"], shuffle=False)
            images, true_labels = next(iter(ldr))
            images = to_device(images)

            initial_net.eval()
            trainer.model.eval()
            with torch.no_grad():
                output_local = initial_net(images).detach()
                output_server = trainer.model(images).detach()
            local_logits = Softmax(output_local / cfg["temp"])
            server_logits = Softmax(output_server / cfg["temp"])
            est_labels, est_idx, est_var, est_ratio = get_label_VAT(
                local_logits, server_logits, cfg["thre"], cfg["comp"])
            total_est_labels += len(est_labels)

            if len(est_labels) != 0:
                ldr_rand = DataLoader(Subset(unsupdataset_rand, indices=data_idx),
                                      batch_size=cfg["bs"], shuffle=False)
                rand_images, _ = next(iter(ldr_rand))
                rand_images = to_device(rand_images)
                est_labels = to_device(est_labels)

                net.train()
                output = net(rand_images[est_idx]) if cfg.get("uda", 0) == 1 \
                    else net(images[est_idx])
                output_norand = net(images[est_idx])

                unsup_loss = loss_func(output, est_labels)
                kl_point = KL_pointLoss(LogSoftmax(output_norand / cfg["temp"]),
                                        LogSoftmax(output_server[est_idx] / cfg["temp"]))
                consist_loss = torch.tensor(0.0, device=output.device)
                consist_tmp = 0
                for i in range(len(est_var)):
                    if torch.argmax(local_logits[est_idx[i]]) == \
                            torch.argmax(server_logits[est_idx[i]]):
                        consist_loss = consist_loss + (kl_point[i] * est_var[i]).sum()
                        consist_tmp += 1
                if consist_tmp:
                    consist_loss = consist_loss / consist_tmp

                reg_loss = torch.tensor(0.0, device=output.device)
                initial_net.eval()
                for p, prev in zip(net.parameters(), initial_net.parameters()):
                    reg_loss = reg_loss + MSELoss(p, prev)

                total = (cfg["unsup_lamb"] * unsup_loss
                         + cfg["vat_consis"] * consist_loss
                         + cfg["l2_lambda"] * reg_loss)
                optimizer.zero_grad()
                total.backward()
                optimizer.step()

    return total_est_labels, sum_train_loss / cfg["ensize"], net.state_dict()

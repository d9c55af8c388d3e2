"""Checkpoint load / resume helpers (reference timm/models/helpers.py:8-93):
`module.` prefix strip, EMA state selection, tolerant resume."""

import logging
import os
from collections import OrderedDict

import torch


def load_state_dict(checkpoint_path, use_ema=False):
    if checkpoint_path and os.path.isfile(checkpoint_path):
        checkpoint = torch.load(checkpoint_path, map_location='cpu',
                                weights_only=False)
        state_dict_key = 'state_dict'
        if isinstance(checkpoint, dict):
            if use_ema and 'state_dict_ema' in checkpoint:
                state_dict_key = 'state_dict_ema'
        if state_dict_key and state_dict_key in checkpoint:
            new_state_dict = OrderedDict()
            for k, v in checkpoint[state_dict_key].items():
                name = k[7:] if k.startswith('module') else k
                new_state_dict[name] = v
            state_dict = new_state_dict
        else:
            state_dict = checkpoint
        logging.info("Loaded %s from checkpoint '%s'", state_dict_key,
                     checkpoint_path)
        return state_dict
    logging.error("No checkpoint found at '%s'", checkpoint_path)
    raise FileNotFoundError()


def load_checkpoint(model, checkpoint_path, use_ema=False, strict=True):
    state_dict = load_state_dict(checkpoint_path, use_ema)
    model.load_state_dict(state_dict, strict=strict)


def resume_checkpoint(model, checkpoint_path):
    other_state = {}
    resume_epoch = None
    if os.path.isfile(checkpoint_path):
        checkpoint = torch.load(checkpoint_path, map_location='cpu',
                                weights_only=False)
        if isinstance(checkpoint, dict) and 'state_dict' in checkpoint:
            new_state_dict = OrderedDict()
            for k, v in checkpoint['state_dict'].items():
                name = k[7:] if k.startswith('module') else k
                new_state_dict[name] = v
            model.load_state_dict(new_state_dict)
            if 'optimizer' in checkpoint:
                other_state['optimizer'] = checkpoint['optimizer']
            if 'amp' in checkpoint:
                other_state['amp'] = checkpoint['amp']
            if 'epoch' in checkpoint:
                resume_epoch = checkpoint['epoch']
                if 'version' in checkpoint and checkpoint['version'] > 1:
                    resume_epoch += 1
            logging.info("Loaded checkpoint '%s' (epoch %s)", checkpoint_path,
                         resume_epoch)
        else:
            model.load_state_dict(checkpoint)
            logging.info("Loaded checkpoint '%s'", checkpoint_path)
        return other_state, resume_epoch
    logging.error("No checkpoint found at '%s'", checkpoint_path)
    raise FileNotFoundError()

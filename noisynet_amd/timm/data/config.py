"""resolve_data_config (reference timm/data/config.py:5-78)."""

import logging

DEFAULT_CROP_PCT = 0.875
IMAGENET_DEFAULT_MEAN = (0., 0., 0.)       # reference normalizes /255 only
IMAGENET_DEFAULT_STD = (255., 255., 255.)


def resolve_data_config(args, default_cfg=None, model=None, verbose=True):
    default_cfg = default_cfg or {}
    if not default_cfg and model is not None and hasattr(model, 'default_cfg'):
        default_cfg = model.default_cfg

    # input size
    in_chans = 3
    if args.get('chk_loc', None):
        pass
    input_size = (in_chans, 224, 224)
    if args.get('img_size') is not None:
        img_size = args['img_size']
        input_size = (in_chans, img_size, img_size)
    elif 'input_size' in default_cfg:
        input_size = default_cfg['input_size']

    data_config = {
        'input_size': input_size,
        'interpolation': args.get('interpolation') or
        default_cfg.get('interpolation', 'bicubic'),
        'mean': args.get('mean') or default_cfg.get('mean', IMAGENET_DEFAULT_MEAN),
        'std': args.get('std') or default_cfg.get('std', IMAGENET_DEFAULT_STD),
        'crop_pct': args.get('crop_pct') or default_cfg.get('crop_pct', DEFAULT_CROP_PCT),
    }
    if verbose:
        logging.info('Data processing configuration: %s', data_config)
    return data_config

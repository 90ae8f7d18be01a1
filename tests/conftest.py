import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: needs a ROCm GPU (run on an MI355X box)')


@pytest.fixture
def base_cfg():
    from heterofl_amd.config import default_config
    cfg = default_config()
    cfg['device'] = 'cpu'
    cfg['engine'] = 'sequential'
    cfg['metric_name'] = {
        'train': {'Local': ['Local-Loss', 'Local-Accuracy']},
        'test': {'Local': ['Local-Loss', 'Local-Accuracy'],
                 'Global': ['Global-Loss', 'Global-Accuracy']}}
    return cfg


def make_cfg(base_cfg, control_name, data_name='MNIST', model_name='conv',
             classes_size=10, num_tokens=64):
    from heterofl_amd.control import process_control, CONTROL_FIELDS
    cfg = dict(base_cfg)
    cfg['control'] = dict(zip(CONTROL_FIELDS, control_name.split('_')))
    cfg['control_name'] = control_name
    cfg['data_name'] = data_name
    cfg['model_name'] = model_name
    process_control(cfg)
    cfg['classes_size'] = classes_size
    if model_name == 'transformer':
        cfg['num_tokens'] = num_tokens
    return cfg

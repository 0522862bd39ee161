import numpy as np
import pytest
import torch

from ding.torch_utils import (
    MLP, fc_block, conv2d_block, one_hot, NoisyLinearLayer, ResBlock, ResFCBlock, LSTM, PytorchLSTM, get_lstm,
    sequence_mask, Transformer, GTrXL, ScatterConnection, PopArt, to_device, to_tensor, to_ndarray, to_list,
    Adam, RMSprop, PCGrad, CheckpointHelper, CountVar, LabelSmoothCELoss, ContrastiveLoss, cov, fold_batch,
    unfold_batch,
)


def test_mlp_and_blocks():
    net = MLP(4, 32, 8, layer_num=3, activation='relu', norm_type='LN')
    x = torch.randn(7, 4)
    assert net(x).shape == (7, 8)
    cb = conv2d_block(3, 16, 3, 1, 1, activation='relu', norm_type='BN')
    assert cb(torch.randn(2, 3, 8, 8)).shape == (2, 16, 8, 8)
    oh = one_hot(torch.tensor([0, 2, -1]), 3)
    assert oh.shape == (3, 3) and oh[2].sum() == 0 and oh[1, 2] == 1


def test_noisy_linear():
    layer = NoisyLinearLayer(8, 4)
    x = torch.randn(5, 8)
    layer.train()
    y1 = layer(x)
    layer.reset_noise()
    y2 = layer(x)
    assert not torch.allclose(y1, y2)
    layer.eval()
    assert torch.allclose(layer(x), layer(x))


def test_res_blocks():
    assert ResBlock(8, norm_type='BN')(torch.randn(2, 8, 6, 6)).shape == (2, 8, 6, 6)
    assert ResBlock(8, norm_type='BN', res_type='downsample')(torch.randn(2, 8, 6, 6)).shape == (2, 8, 3, 3)
    assert ResFCBlock(16)(torch.randn(3, 16)).shape == (3, 16)


@pytest.mark.parametrize('lstm_type', ['normal', 'pytorch', 'gru'])
def test_lstm_variants(lstm_type):
    T, B, C, H = 4, 3, 8, 16
    net = get_lstm(lstm_type, C, H, num_layers=2)
    x = torch.randn(T, B, C)
    out, state = net(x, None, list_next_state=True)
    assert out.shape == (T, B, H)
    assert len(state) == B and 'h' in state[0]
    out2, state2 = net(x, state, list_next_state=False)
    assert out2.shape == (T, B, H)
    # grad flows
    out2.mean().backward()


def test_lstm_statefulness():
    net = LSTM(4, 8, 1)
    x = torch.randn(2, 1, 4)
    _, s1 = net(x, None, list_next_state=False)
    out_a, _ = net(x, s1, list_next_state=False)
    out_b, _ = net(x, None, list_next_state=False)
    assert not torch.allclose(out_a, out_b)


def test_sequence_mask():
    m = sequence_mask(torch.tensor([1, 3]), max_len=4)
    assert m.tolist() == [[True, False, False, False], [True, True, True, False]]


def test_transformer():
    net = Transformer(input_dim=16, head_dim=8, hidden_dim=32, output_dim=32, head_num=2, mlp_num=2, layer_num=2)
    x = torch.randn(2, 5, 16)
    mask = torch.ones(2, 5, dtype=torch.bool)
    assert net(x, mask).shape == (2, 5, 32)


def test_gtrxl_memory():
    net = GTrXL(input_dim=12, head_dim=8, embedding_dim=16, head_num=2, mlp_num=2, layer_num=2, memory_len=6)
    x = torch.randn(4, 3, 12)
    out = net(x)
    assert out['logit'].shape == (4, 3, 16)
    assert out['memory'].shape == (3, 6, 3, 16)  # [L+1, mem, B, C]
    out2 = net(torch.randn(4, 3, 12))
    assert not torch.allclose(out['memory'], out2['memory'])


def test_scatter_connection():
    sc = ScatterConnection('add')
    B, M, N, H, W = 2, 4, 3, 5, 6
    x = torch.randn(B, M, N)
    loc = torch.stack([torch.randint(0, H, (B, M)), torch.randint(0, W, (B, M))], dim=-1)
    out = sc(x, (H, W), loc)
    assert out.shape == (B, N, H, W)
    assert torch.allclose(out.sum(), x.sum(), atol=1e-5)
    sc2 = ScatterConnection('cover')
    assert sc2(x, (H, W), loc).shape == (B, N, H, W)


def test_popart():
    head = PopArt(8, 1)
    x = torch.randn(16, 8)
    out = head(x)
    before = head(x)['unnormalized_pred']
    head.update_parameters(torch.randn(16, 1) * 10 + 5)
    after = head(x)['unnormalized_pred']
    assert torch.allclose(before, after, atol=1e-3)  # output-preserving


def test_data_helpers():
    d = {'a': np.ones((2, 2)), 'b': [1, 2], 'c': 1.5}
    t = to_tensor(d)
    assert isinstance(t['a'], torch.Tensor) and t['a'].dtype == torch.float32
    n = to_ndarray(t)
    assert isinstance(n['a'], np.ndarray)
    l = to_list(t)
    assert l['b'] == [1.0, 2.0]
    assert to_device(t, 'cpu')['a'].device.type == 'cpu'


def test_optimizers():
    net = torch.nn.Linear(4, 2)
    opt = Adam(net.parameters(), lr=1e-3, grad_clip_type='clip_norm', clip_value=0.5)
    loss = net(torch.randn(8, 4)).pow(2).mean() * 1000
    loss.backward()
    opt.step()
    from ding.torch_utils import calculate_grad_norm
    net2 = torch.nn.Linear(4, 2)
    opt2 = RMSprop(net2.parameters(), lr=1e-3, grad_clip_type='ignore_norm', ignore_value=1e-9)
    loss2 = net2(torch.randn(8, 4)).pow(2).mean() * 1000
    loss2.backward()
    w_before = net2.weight.clone()
    opt2.step()
    assert torch.allclose(net2.weight, w_before)  # grads zeroed by ignore


def test_pcgrad():
    net = torch.nn.Linear(4, 2)
    opt = PCGrad(torch.optim.SGD(net.parameters(), lr=0.1))
    x = torch.randn(8, 4)
    losses = [net(x)[:, 0].mean(), -net(x)[:, 0].mean() + net(x)[:, 1].mean()]
    opt.pc_backward(losses)
    opt.step()


def test_checkpoint_helper(tmp_path):
    net = torch.nn.Linear(3, 3)
    helper = CheckpointHelper()
    it = CountVar(7)
    p = str(tmp_path / "ckpt.pth.tar")
    helper.save(p, net, optimizer=torch.optim.Adam(net.parameters()), last_iter=it)
    net2 = torch.nn.Linear(3, 3)
    it2 = CountVar(0)
    helper.load(p, net2, last_iter=it2)
    assert it2.val == 7
    assert torch.equal(net.weight, net2.weight)


def test_losses_and_math():
    ce = LabelSmoothCELoss(0.1)
    loss = ce(torch.randn(4, 5), torch.tensor([0, 1, 2, 3]))
    assert loss.shape == ()
    cl = ContrastiveLoss(8, 8)
    assert cl(torch.randn(6, 8), torch.randn(6, 8)).shape == ()
    x = torch.randn(10, 3)
    c = cov(x)
    assert np.allclose(c.numpy(), np.cov(x.numpy(), rowvar=False), atol=1e-5)
    folded, dims = fold_batch(torch.randn(4, 3, 7))
    assert folded.shape == (12, 7)
    assert unfold_batch(folded, dims).shape == (4, 3, 7)

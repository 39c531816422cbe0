"""Model zoo forward-shape tests (CPU)."""
import pytest
import torch

from mgwfbp_amd import models

CIFAR = ['resnet20', 'resnet56', 'resnet110', 'resnet_mod20',
         'preresnet20', 'preresnet110', 'resnext29_8_64',
         'densenet100_12', 'caffe_cifar', 'vgg16', 'vgg19']
IMAGENET = ['resnet18', 'resnet50', 'alexnet', 'googlenet', 'vgg16i']


@pytest.mark.parametrize('name', CIFAR)
def test_cifar_models(name):
    net = getattr(models, name)()
    y = net(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)


@pytest.mark.parametrize('name', IMAGENET)
def test_imagenet_models(name):
    net = getattr(models, name)()
    y = net(torch.randn(1, 3, 224, 224))
    assert y.shape == (1, 1000)


def test_resnet50_param_count():
    # canonical ResNet-50 = 25,557,032 parameters
    net = models.resnet50()
    assert sum(p.numel() for p in net.parameters()) == 25557032


def test_inception_models():
    assert models.inceptionv3()(torch.randn(1, 3, 299, 299)).shape == \
        (1, 1000)
    assert models.inceptionv4()(torch.randn(1, 3, 299, 299)).shape == \
        (1, 1000)


def test_small_models():
    assert models.LeNet()(torch.randn(2, 3, 32, 32)).shape == (2, 10)
    assert models.MnistNet()(torch.randn(2, 1, 28, 28)).shape == (2, 10)
    assert models.FCN5Net()(torch.randn(2, 1, 28, 28)).shape == (2, 10)
    assert models.LinearRegression()(torch.randn(2, 1, 28, 28)).shape == (2, 10)


def test_ptb_lstm():
    m = models.lstm(vocab_size=1000, batch_size=2, embedding_dim=64,
                    hidden_dim=64)
    h = m.init_hidden()
    out, h2 = m(torch.randint(0, 1000, (2, 35)), h)
    assert out.shape == (2, 35, 1000)
    h3 = models.repackage_hidden(h2)
    assert not h3[0].requires_grad


def test_deepspeech_lengths():
    net, ext = models.LSTMAN4(rnn_hidden_size=64, nb_layers=2)
    x = torch.randn(2, 1, 161, 120)
    lens = torch.tensor([120, 80])
    out, olens = net(x, lens)
    assert out.shape[1] == 2 and out.shape[2] == 29
    assert olens[0] == out.shape[0]          # T' of the longest
    assert olens[1] < olens[0]


def test_greedy_decoder_collapses_repeats_and_blanks():
    dec = models.GreedyDecoder()
    # logits favoring _ A A _ B  -> "AB"
    T, C = 5, 29
    logits = torch.full((1, T, C), -10.0)
    # LABELS = "_'ABC..." — blank=0, repeats collapse: _ ' ' _ A -> "'A"
    seq = [0, 1, 1, 0, 2]
    for t, c in enumerate(seq):
        logits[0, t, c] = 10.0
    out = dec.decode(logits)
    assert out[0] == "'A"


def test_wer():
    dec = models.GreedyDecoder()
    assert dec.wer('a b c', 'a b c') == 0
    assert dec.wer('a x c', 'a b c') == pytest.approx(1 / 3)

"""The examples train (parity with the reference's examples/mnist, iris)."""
import runpy
import sys


def test_iris_example():
    sys.path.insert(0, "examples")
    from examples.iris import main
    assert main(steps=120) > 0.9


def test_mnist_example():
    from examples.mnist_ffnn import main
    acc = main(epochs=2)
    assert acc > 0.3   # well above 10% chance on 10 classes

"""The reference's small e2e workload families (test/mnist/*.yaml,
test/tensorflow/t1.yaml LSTM, test/cifar10/): a compact CNN and an
LSTM classifier. They exist for workload parity — fractional pods in
the reference's own tests are mostly these small jobs, which exercise
the bursty/idle-release side of the token scheduler rather than the
saturating side ResNet50 covers."""
import torch
import torch.nn as nn


class MnistCNN(nn.Module):
    """Conv-Conv-FC digit classifier (28x28x1 inputs)."""

    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(1, 32, 3, padding=1), nn.ReLU(inplace=True),
            nn.MaxPool2d(2),
            nn.Conv2d(32, 64, 3, padding=1), nn.ReLU(inplace=True),
            nn.MaxPool2d(2),
        )
        self.classifier = nn.Sequential(
            nn.Flatten(), nn.Linear(64 * 7 * 7, 128),
            nn.ReLU(inplace=True), nn.Linear(128, num_classes),
        )

    def forward(self, x):
        return self.classifier(self.features(x))


class LSTMClassifier(nn.Module):
    """Sequence classifier (the reference's t1.yaml LSTM workload
    shape: batch of sequences -> hidden state -> logits)."""

    def __init__(self, vocab: int = 1000, embed: int = 128,
                 hidden: int = 256, num_classes: int = 10,
                 layers: int = 2):
        super().__init__()
        self.embedding = nn.Embedding(vocab, embed)
        self.lstm = nn.LSTM(embed, hidden, num_layers=layers,
                            batch_first=True)
        self.head = nn.Linear(hidden, num_classes)

    def forward(self, tokens):
        x = self.embedding(tokens)
        out, _ = self.lstm(x)
        return self.head(out[:, -1])


def mnist_cnn(num_classes: int = 10):
    return MnistCNN(num_classes)


def lstm(num_classes: int = 10):
    return LSTMClassifier(num_classes=num_classes)


def synthetic_batch(model: nn.Module, batch: int = 32,
                    device: str = "cpu"):
    """Matching (input, target) synthetic batch for either family."""
    if isinstance(model, LSTMClassifier):
        x = torch.randint(0, model.embedding.num_embeddings,
                          (batch, 64), device=device)
        y = torch.randint(0, model.head.out_features, (batch,),
                          device=device)
    else:
        x = torch.randn(batch, 1, 28, 28, device=device)
        y = torch.randint(0, 10, (batch,), device=device)
    return x, y

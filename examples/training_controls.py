"""Training controls: EarlyStopping callbacks, validation_split and the
engine's val_loss history — the keras-config surface on the MI355X
engine.

Run: python examples/training_controls.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from gordo_amd.machine.model.models import KerasAutoEncoder


def main():
    rng = np.random.default_rng(0)
    X = rng.random((512, 16))

    model = KerasAutoEncoder(
        kind="feedforward_hourglass",
        epochs=100,                  # ceiling; EarlyStopping decides
        batch_size=64,
        validation_split=0.2,        # keras semantics: last 20% held out
        callbacks=[{
            "tensorflow.keras.callbacks.EarlyStopping": {
                "monitor": "val_loss",
                "patience": 3,
                "min_delta": 1e-3,
            }
        }],
    )
    model.fit(X)
    hist = model.get_metadata()["history"]
    print(f"stopped after {len(hist['loss'])} epochs (ceiling was 100)")
    print(f"final loss {hist['loss'][-1]:.5f}  "
          f"val_loss {hist['val_loss'][-1]:.5f}")
    assert len(hist["loss"]) < 100
    assert np.isfinite(hist["val_loss"]).all()


if __name__ == "__main__":
    main()

"""Importable serving app for the MNIST-shape MLP (generalized kernels).

`unionml-amd serve unionml_amd.models.mnist_serve:app --model-path M` —
usable with `--workers N` for multi-process serving (each worker loads
the artifact from $UNIONML_MODEL_PATH and, on an MI355X, captures its
own bucketed inference hipGraphs over the generalized predict kernel
at startup).
"""

import torch
from fastapi import FastAPI

from unionml_amd.models.mnist import model

app = FastAPI()
model.serve(app, batch=torch.cuda.is_available())

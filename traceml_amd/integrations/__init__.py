"""Framework integrations: HuggingFace Trainer, PyTorch Lightning, Ray
Train, Accelerate. Each adapter declares the step-time streams it owes
(see _capability) and is covered by the stream-conformance test gate."""

"""Reference-layout alias package (Theano-MPI wrapped a Keras Wide-ResNet
under models/keras_model_zoo/ to prove the model contract is
framework-agnostic — SURVEY.md §2.4)."""

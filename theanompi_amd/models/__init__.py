"""Model zoo + the model contract.

The model duck-type contract (Theano-MPI README / SURVEY.md §1 L4->L2 —
"the single most important API to preserve"): a model class takes a
``config`` dict (rank/size/device/verbose ...) and exposes

  attributes : params (list of trainable tensors), data (data object),
               batch_size, n_epochs, current_info fields
  methods    : compile_iter_fns(sync_type), train_iter(count, recorder),
               val_iter(count, recorder), adjust_hyperp(epoch),
               load(path, epoch), save(path, epoch), reset_iter(mode)

Any object satisfying it trains under any rule (BSP/EASGD/GOSGD) — the
reference proved the point by wrapping Keras and Lasagne models
(SURVEY.md §2.4); here the zoo models share a common base class
(models/base_model.py) but the workers only touch the contract.
"""

"""Reference-layout alias package (Theano-MPI kept Lasagne-recipe models
under models/lasagne_model_zoo/ — SURVEY.md §2.4).  Here the same
architectures are native citizens of the unified layer library; these
modules re-export them so reference session scripts that import
theanompi.models.lasagne_model_zoo.* keep working."""

// Runtime bindings: planner / scheduler / executor / MPI / state / snapshot.
// Filled in as each subsystem lands.
#include <pybind11/pybind11.h>

namespace py = pybind11;

void initRuntimeBindings(py::module_& m)
{
    (void)m;
}

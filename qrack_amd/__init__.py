"""qrack_amd — MI355X-native quantum computer simulator.

A from-scratch framework with the capabilities of unitaryfoundation/qrack
(QInterface API, layer stack, save/load formats), built for AMD MI355X
(gfx950): hand-written HIP/CDNA4 kernels for the state-vector hot path and
torch.distributed (RCCL over xGMI) for multi-GPU paging.

Reference parity map (see SURVEY.md at repo root): the Python-visible
simulator class corresponds to the reference's pinvoke simulator handle
(/root/reference/src/pinvoke_api.cpp), and layer names to the
QInterfaceEngine enum (/root/reference/include/qinterface.hpp:37-132).
"""

from qrack_amd._qrack import (  # noqa: F401
    QInterfaceF,
    QInterfaceD,
    QCircuitF,
    QCircuitD,
    QNeuronF,
    QNeuronD,
    create,
    create_d,
    hip_device_count,
    profile_report,
    profile_reset,
    save_stabilizer_F,
    save_stabilizer_D,
    load_stabilizer_F,
    load_stabilizer_D,
    lossy_save_F,
    lossy_save_D,
    lossy_load_F,
    lossy_load_D,
    __version__,
)

Pauli_I, Pauli_X, Pauli_Z, Pauli_Y = 0, 1, 2, 3


def create_simulator(
    qubits,
    precision="fp32",
    layers=None,
    engine="auto",
    init_perm=0,
    seed=-1,
    device_id=-1,
    pages_per_device=1,
    devices=None,
):
    """Create a simulator stack.

    engine="auto" picks "hip" when a GPU is visible, else "cpu".
    layers, when given, is the explicit outer-to-inner layer list
    (e.g. ["pager", "hip"]).
    """
    if layers == "optimal":
        # the canonical arranged stack (reference: CreateArrangedLayersFull,
        # qfactory.hpp:266-314): Schmidt decomposition over tableau-hybrid
        # over the width-switched CPU/GPU engine
        layers = ["qunit", "stabilizer_hybrid", "hybrid"]
    if layers is None:
        if engine == "auto":
            engine = "hip" if hip_device_count() > 0 else "cpu"
        layers = [engine]
    fn = create if precision == "fp32" else create_d
    return fn(
        qubits,
        layers=list(layers),
        init_perm=init_perm,
        seed=seed,
        device_id=device_id,
        pages_per_device=pages_per_device,
        devices=list(devices) if devices else [],
    )

from .sh import (
    get_spherical_harmonics,
    get_spherical_harmonics_element,
    clear_spherical_harmonics_cache,
    sh_packed_from_angles,
    sh_packed_from_cartesian,
)
from .wigner import rot, rot_y, rot_z, compose, irr_repr, spherical_harmonics, x_to_alpha_beta, wigner_d
from .basis import basis_transformation_Q_J, get_basis, get_basis_packed, get_R_tensor

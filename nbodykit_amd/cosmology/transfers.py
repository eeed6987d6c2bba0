"""
Eisenstein & Hu (1998) matter transfer functions (fitting formulas from
the published paper, "Baryonic Features in the Matter Transfer Function",
ApJ 496, 605).  Restates ``nbodykit/cosmology/power/transfers.py:73-255``:
normalized to unity as k -> 0 at z = 0, with redshift scaling through
``Cosmology.scale_independent_growth_factor``.
"""
import numpy


class EisensteinHu(object):
    """EH98 transfer with BAO wiggles (paper eqs. 2-24; reference
    transfers.py:73-182)."""

    def __init__(self, cosmo, redshift):
        self.cosmo = cosmo
        self.redshift = redshift

        Obh2 = cosmo.Omega0_b * cosmo.h ** 2
        Omh2 = cosmo.Omega0_m * cosmo.h ** 2
        fb = cosmo.Omega0_b / cosmo.Omega0_m
        theta = cosmo.Tcmb0 / 2.7
        self.Obh2, self.Omh2, self.f_baryon, self.theta_cmb = Obh2, Omh2, fb, theta

        # equality epoch (EH98 eqs. 2-3); z_eq here is 1+z_eq
        self.z_eq = 2.5e4 * Omh2 * theta ** -4
        self.k_eq = 0.0746 * Omh2 * theta ** -2          # 1/Mpc

        # drag epoch (eq. 4)
        b1 = 0.313 * Omh2 ** -0.419 * (1 + 0.607 * Omh2 ** 0.674)
        b2 = 0.238 * Omh2 ** 0.223
        self.z_drag = (1291 * Omh2 ** 0.251 / (1. + 0.659 * Omh2 ** 0.828)
                       * (1. + b1 * Obh2 ** b2))

        # baryon-to-photon momentum ratio (eq. 5) and sound horizon (eq. 6)
        self.r_drag = 31.5 * Obh2 * theta ** -4 * (1000. / (1 + self.z_drag))
        self.r_eq = 31.5 * Obh2 * theta ** -4 * (1000. / self.z_eq)
        self.sound_horizon = (2. / (3. * self.k_eq)
                              * numpy.sqrt(6. / self.r_eq)
                              * numpy.log((numpy.sqrt(1 + self.r_drag)
                                           + numpy.sqrt(self.r_drag + self.r_eq))
                                          / (1 + numpy.sqrt(self.r_eq))))

        # Silk damping scale (eq. 7)
        self.k_silk = (1.6 * Obh2 ** 0.52 * Omh2 ** 0.73
                       * (1 + (10.4 * Omh2) ** -0.95))

        # CDM suppression alpha_c (eq. 11)
        a1 = (46.9 * Omh2) ** 0.670 * (1 + (32.1 * Omh2) ** -0.532)
        a2 = (12.0 * Omh2) ** 0.424 * (1 + (45.0 * Omh2) ** -0.582)
        self.alpha_c = a1 ** -fb * a2 ** (-fb ** 3)

        # CDM shift beta_c (eq. 12)
        bc1 = 0.944 / (1 + (458 * Omh2) ** -0.708)
        bc2 = 0.395 * Omh2 ** -0.0266
        self.beta_c = 1. / (1 + bc1 * ((1 - fb) ** bc2) - 1)

        # baryon envelope alpha_b (eqs. 14-15)
        y = self.z_eq / (1 + self.z_drag)
        G = y * (-6. * numpy.sqrt(1 + y)
                 + (2. + 3. * y) * numpy.log((numpy.sqrt(1 + y) + 1)
                                             / (numpy.sqrt(1 + y) - 1)))
        self.alpha_b = (2.07 * self.k_eq * self.sound_horizon
                        * (1 + self.r_drag) ** -0.75 * G)

        # baryon oscillation node shift (eqs. 23-24)
        self.beta_node = 8.41 * Omh2 ** 0.435
        self.beta_b = (0.5 + fb
                       + (3. - 2. * fb) * numpy.sqrt((17.2 * Omh2) ** 2 + 1))

    def __call__(self, k):
        """T(k) for k in h/Mpc, normalized to 1 at k->0, scaled by D1(z)."""
        if numpy.isscalar(k) and k == 0.:
            return 1.0

        k = numpy.asarray(k)
        valid = k > 0.

        kMpc = k[valid] * self.cosmo.h     # 1/Mpc
        q = kMpc / (13.41 * self.k_eq)
        ks = kMpc * self.sound_horizon

        # CDM piece (eqs. 17-20)
        ln_beta = numpy.log(numpy.e + 1.8 * self.beta_c * q)
        ln_nobeta = numpy.log(numpy.e + 1.8 * q)
        C_alpha = 14.2 / self.alpha_c + 386. / (1 + 69.9 * q ** 1.08)
        C_noalpha = 14.2 + 386. / (1 + 69.9 * q ** 1.08)

        def T0(a, b):
            return a / (a + b * q ** 2)

        f = 1. / (1. + (ks / 5.4) ** 4)
        T_c = f * T0(ln_beta, C_noalpha) + (1 - f) * T0(ln_beta, C_alpha)

        # baryon piece (eqs. 21-24)
        s_tilde = self.sound_horizon * (1 + (self.beta_node / ks) ** 3) ** (-1. / 3.)
        T_b = (T0(ln_nobeta, C_noalpha) / (1 + (ks / 5.2) ** 2)
               + self.alpha_b / (1 + (self.beta_b / ks) ** 3)
               * numpy.exp(-(kMpc / self.k_silk) ** 1.4))
        T_b = numpy.sinc(kMpc * s_tilde / numpy.pi) * T_b

        T = numpy.ones(valid.shape)
        T[valid] = self.f_baryon * T_b + (1 - self.f_baryon) * T_c
        return T * self.cosmo.scale_independent_growth_factor(self.redshift)


class NoWiggleEisensteinHu(object):
    """EH98 zero-baryon ("no-wiggle") shape fit (paper eqs. 28-31;
    reference transfers.py:184-255)."""

    def __init__(self, cosmo, redshift):
        self.cosmo = cosmo
        self.redshift = redshift

        Obh2 = cosmo.Omega0_b * cosmo.h ** 2
        Omh2 = cosmo.Omega0_m * cosmo.h ** 2
        fb = cosmo.Omega0_b / cosmo.Omega0_m
        theta = cosmo.Tcmb0 / 2.7
        self.Obh2, self.Omh2, self.f_baryon, self.theta_cmb = Obh2, Omh2, fb, theta

        self.k_eq = 0.0746 * Omh2 * theta ** -2          # 1/Mpc
        # approximate sound horizon (eq. 26), in Mpc/h
        self.sound_horizon = (cosmo.h * 44.5 * numpy.log(9.83 / Omh2)
                              / numpy.sqrt(1 + 10 * Obh2 ** 0.75))
        # shape-parameter suppression (eq. 31)
        self.alpha_gamma = (1 - 0.328 * numpy.log(431 * Omh2) * fb
                            + 0.38 * numpy.log(22.3 * Omh2) * fb ** 2)

    def __call__(self, k):
        if numpy.isscalar(k) and k == 0.:
            return 1.0

        k = numpy.asarray(k)
        valid = k > 0.

        kMpc = k[valid] * self.cosmo.h      # 1/Mpc
        ks = kMpc * self.sound_horizon / self.cosmo.h
        q = kMpc / (13.41 * self.k_eq)

        gamma_eff = self.Omh2 * (self.alpha_gamma
                                 + (1 - self.alpha_gamma) / (1 + (0.43 * ks) ** 4))
        q_eff = q * self.Omh2 / gamma_eff
        L0 = numpy.log(2 * numpy.e + 1.8 * q_eff)
        C0 = 14.2 + 731.0 / (1 + 62.5 * q_eff)

        T = numpy.ones(valid.shape)
        T[valid] = L0 / (L0 + C0 * q_eff ** 2)
        return T * self.cosmo.scale_independent_growth_factor(self.redshift)

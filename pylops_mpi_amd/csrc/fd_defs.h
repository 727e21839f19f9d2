// fd_defs.h — shared finite-difference stencil tables, halo accessor and
// edge fixups used by both the distributed (pam.hip) and serial
// (fdserial.hip) kernels.  See pam.hip header comment for design notes.
#ifndef PAM_FD_DEFS_H
#define PAM_FD_DEFS_H

#include <hip/hip_runtime.h>
#include <stdint.h>

template <typename T> struct VecW;                // 16-B vector width per T
template <> struct VecW<double> { static constexpr int value = 2; };
template <> struct VecW<float> { static constexpr int value = 4; };

template <typename T, int V> struct VecT;
template <> struct VecT<double, 2> { using type = double2; };
template <> struct VecT<double, 4> { using type = double4; };
template <> struct VecT<float, 2> { using type = float2; };
template <> struct VecT<float, 4> { using type = float4; };

template <typename T, int V>
__device__ __forceinline__ void loadv(const T* __restrict__ p, T* v) {
  if constexpr (V == 1) {
    v[0] = *p;
  } else if constexpr (sizeof(T) == 8 && V == 4) {
    // two 16-B loads (32 B/lane; no native 32-B vector load)
    double2 a = *reinterpret_cast<const double2*>(p);
    double2 b = *reinterpret_cast<const double2*>(p + 2);
    v[0] = a.x;
    v[1] = a.y;
    v[2] = b.x;
    v[3] = b.y;
  } else {
    using VT = typename VecT<T, V>::type;
    VT t = *reinterpret_cast<const VT*>(p);
    v[0] = t.x;
    v[1] = t.y;
    if constexpr (V == 4) {
      v[2] = t.z;
      v[3] = t.w;
    }
  }
}

// nontemporal-load variant (TH_NT): the rolling stencil reads every
// input element exactly once, so L2 allocation on its loads is pure
// overhead — A/B'd via PAM_FD_NTL (r02).
template <typename T, int V>
__device__ __forceinline__ void loadv_nt(const T* __restrict__ p, T* v) {
  if constexpr (V == 1) {
    v[0] = __builtin_nontemporal_load(p);
  } else if constexpr (sizeof(T) == 8 && V == 4) {
    v[0] = __builtin_nontemporal_load(p);
    v[1] = __builtin_nontemporal_load(p + 1);
    v[2] = __builtin_nontemporal_load(p + 2);
    v[3] = __builtin_nontemporal_load(p + 3);
  } else {
#pragma unroll
    for (int k = 0; k < V; ++k) v[k] = __builtin_nontemporal_load(p + k);
  }
}

template <typename T, int V>
__device__ __forceinline__ void storev(T* __restrict__ p, const T* v) {
  if constexpr (V == 1) {
    *p = v[0];
  } else if constexpr (sizeof(T) == 8 && V == 4) {
    double2 a, b;
    a.x = v[0];
    a.y = v[1];
    b.x = v[2];
    b.y = v[3];
    *reinterpret_cast<double2*>(p) = a;
    *reinterpret_cast<double2*>(p + 2) = b;
  } else {
    using VT = typename VecT<T, V>::type;
    VT t;
    t.x = v[0];
    t.y = v[1];
    if constexpr (V == 4) {
      t.z = v[2];
      t.w = v[3];
    }
    *reinterpret_cast<VT*>(p) = t;
  }
}

// non-temporal (nt cache hint) store for the streaming stencil output.
// Measured (r01): +9% in the standalone probe (scripts/probe_nt_store.hip,
// weak launch geometry) but NO change on the production kernel, which is
// already at the read+write mix ceiling — kept behind PAM_FD_NT=1 as a
// documented negative; nt LOADS are -25% (row re-use lives in L2).
typedef double __attribute__((ext_vector_type(2))) ntv_d2;
typedef float __attribute__((ext_vector_type(4))) ntv_f4;
template <typename T> struct NtVec;
template <> struct NtVec<double> { using type = ntv_d2; static constexpr int W = 2; };
template <> struct NtVec<float> { using type = ntv_f4; static constexpr int W = 4; };

template <typename T, int V>
__device__ __forceinline__ void storev_nt(T* __restrict__ p, const T* v) {
  if constexpr (V < NtVec<T>::W) {  // incl. V==1 and PAM_FD_VEC=2 float
#pragma unroll
    for (int k = 0; k < V; ++k) __builtin_nontemporal_store(v[k], p + k);
  } else {
    // 16-B native-vector nt stores (clang ext_vector: the builtin rejects
    // HIP_vector_type)
    using VT = typename NtVec<T>::type;
    constexpr int W = NtVec<T>::W;
#pragma unroll
    for (int c = 0; c < V / W; ++c) {
      VT t;
#pragma unroll
      for (int k = 0; k < W; ++k) t[k] = v[c * W + k];
      __builtin_nontemporal_store(t, reinterpret_cast<VT*>(p) + c);
    }
  }
}

template <typename T, int V, bool NT>
__device__ __forceinline__ void storev_p(T* __restrict__ p, const T* v) {
  if constexpr (NT)
    storev_nt<T, V>(p, v);
  else
    storev<T, V>(p, v);
}


struct Term {
  int off;       // row offset of the input sample
  double coeff;  // stencil coefficient
  int lo;        // active iff lo <= g <= N-1-hi
  int hi;
};

template <int OP> struct FDDef;
// fd1 forward matvec: y_g = (x_{g+1} - x_g),  g <= N-2      (ref :141-150)
template <> struct FDDef<0> {
  static constexpr int NT = 2, W = 1;
  static constexpr Term TERMS[NT] = {{1, 1.0, 0, 1}, {0, -1.0, 0, 1}};
};
// fd1 forward rmatvec: y_g = x_{g-1}[g>=1] - x_g[g<=N-2]    (ref :153-168)
template <> struct FDDef<1> {
  static constexpr int NT = 2, W = 1;
  static constexpr Term TERMS[NT] = {{0, -1.0, 0, 1}, {-1, 1.0, 1, 0}};
};
// fd1 backward matvec: y_g = (x_g - x_{g-1}), g >= 1        (ref :171-180)
template <> struct FDDef<2> {
  static constexpr int NT = 2, W = 1;
  static constexpr Term TERMS[NT] = {{0, 1.0, 1, 0}, {-1, -1.0, 1, 0}};
};
// fd1 backward rmatvec: y_g = -x_{g+1}[g<=N-2] + x_g[g>=1]  (ref :183-198)
template <> struct FDDef<3> {
  static constexpr int NT = 2, W = 1;
  static constexpr Term TERMS[NT] = {{1, -1.0, 0, 1}, {0, 1.0, 1, 0}};
};
// fd1 centered3 matvec: y_g = 0.5(x_{g+1}-x_{g-1}), 1<=g<=N-2 (ref :201-218)
template <> struct FDDef<4> {
  static constexpr int NT = 2, W = 1;
  static constexpr Term TERMS[NT] = {{1, 0.5, 1, 1}, {-1, -0.5, 1, 1}};
};
// fd1 centered3 rmatvec: y_g = -0.5 x_{g+1}[g<=N-3] + 0.5 x_{g-1}[g>=2]
//                                                            (ref :221-246)
template <> struct FDDef<5> {
  static constexpr int NT = 2, W = 1;
  static constexpr Term TERMS[NT] = {{1, -0.5, 0, 2}, {-1, 0.5, 2, 0}};
};
// fd1 centered5 matvec, 2<=g<=N-3                            (ref :249-273)
template <> struct FDDef<6> {
  static constexpr int NT = 4, W = 2;
  static constexpr Term TERMS[NT] = {{-2, 1.0 / 12, 2, 2},
                                     {-1, -2.0 / 3, 2, 2},
                                     {1, 2.0 / 3, 2, 2},
                                     {2, -1.0 / 12, 2, 2}};
};
// fd1 centered5 rmatvec                                      (ref :276-318)
template <> struct FDDef<7> {
  static constexpr int NT = 4, W = 2;
  static constexpr Term TERMS[NT] = {{2, 1.0 / 12, 0, 4},
                                     {1, -2.0 / 3, 1, 3},
                                     {-1, 2.0 / 3, 3, 1},
                                     {-2, -1.0 / 12, 4, 0}};
};
// fd2 forward matvec: y_g = x_{g+2}-2x_{g+1}+x_g, g<=N-3     (ref fd2 :124-133)
template <> struct FDDef<8> {
  static constexpr int NT = 3, W = 2;
  static constexpr Term TERMS[NT] = {{2, 1.0, 0, 2},
                                     {1, -2.0, 0, 2},
                                     {0, 1.0, 0, 2}};
};
// fd2 forward rmatvec                                        (ref fd2 :135-160)
template <> struct FDDef<9> {
  static constexpr int NT = 3, W = 2;
  static constexpr Term TERMS[NT] = {{0, 1.0, 0, 2},
                                     {-1, -2.0, 1, 1},
                                     {-2, 1.0, 2, 0}};
};
// fd2 backward matvec: y_g = x_g-2x_{g-1}+x_{g-2}, g>=2      (ref fd2 :162-172)
template <> struct FDDef<10> {
  static constexpr int NT = 3, W = 2;
  static constexpr Term TERMS[NT] = {{0, 1.0, 2, 0},
                                     {-1, -2.0, 2, 0},
                                     {-2, 1.0, 2, 0}};
};
// fd2 backward rmatvec                                       (ref fd2 :174-199)
template <> struct FDDef<11> {
  static constexpr int NT = 3, W = 2;
  static constexpr Term TERMS[NT] = {{2, 1.0, 0, 2},
                                     {1, -2.0, 1, 1},
                                     {0, 1.0, 2, 0}};
};
// fd2 centered matvec: y_g = x_{g+1}-2x_g+x_{g-1}, 1<=g<=N-2 (ref fd2 :201-219)
// W=2: the edge branch reads offsets +-2 (ref fd2 :213-217).
template <> struct FDDef<12> {
  static constexpr int NT = 3, W = 2;
  static constexpr Term TERMS[NT] = {{1, 1.0, 1, 1},
                                     {0, -2.0, 1, 1},
                                     {-1, 1.0, 1, 1}};
};
// fd2 centered rmatvec                                       (ref fd2 :221-256)
template <> struct FDDef<13> {
  static constexpr int NT = 3, W = 2;
  static constexpr Term TERMS[NT] = {{1, 1.0, 0, 2},
                                     {0, -2.0, 1, 1},
                                     {-1, 1.0, 2, 0}};
};


template <typename T>
struct Rows {
  const T* __restrict__ x;
  const T* __restrict__ gf;  // [w, m] trailing planes of rank-1 (or null)
  const T* __restrict__ gb;  // [w, m] leading planes of rank+1 (or null)
  int64_t nloc, m;
  int w;
  __device__ __forceinline__ const T* row(int64_t i) const {
    if (i < 0) return gf + (i + w) * m;
    if (i >= nloc) return gb + (i - nloc) * m;
    return x + i * m;
  }
};

// edge fixups (ref FirstDerivative.py:212-216,238-244,265-271,308-316;
// SecondDerivative.py:213-217,246-254).  Offsets are relative to row i and
// stay within the op's halo width.
template <typename T, int OP, int V>
__device__ __forceinline__ void fd_edge(const Rows<T>& R, int64_t i, int64_t j,
                                        int64_t g, int64_t N, T* acc) {
  T u[V], v[V], w_[V];
  if constexpr (OP == 4) {  // c3 matvec: overwrite boundary rows
    if (g == 0) {
      loadv<T, V>(R.row(i + 1) + j, u);
      loadv<T, V>(R.row(i) + j, v);
      for (int k = 0; k < V; ++k) acc[k] = u[k] - v[k];
    } else if (g == N - 1) {
      loadv<T, V>(R.row(i) + j, u);
      loadv<T, V>(R.row(i - 1) + j, v);
      for (int k = 0; k < V; ++k) acc[k] = u[k] - v[k];
    }
  } else if constexpr (OP == 5) {  // c3 rmatvec: additive
    if (g == 0) {
      loadv<T, V>(R.row(i) + j, u);
      for (int k = 0; k < V; ++k) acc[k] -= u[k];
    } else if (g == 1) {
      loadv<T, V>(R.row(i - 1) + j, u);
      for (int k = 0; k < V; ++k) acc[k] += u[k];
    }
    if (g == N - 2) {
      loadv<T, V>(R.row(i + 1) + j, u);
      for (int k = 0; k < V; ++k) acc[k] -= u[k];
    } else if (g == N - 1) {
      loadv<T, V>(R.row(i) + j, u);
      for (int k = 0; k < V; ++k) acc[k] += u[k];
    }
  } else if constexpr (OP == 6) {  // c5 matvec: overwrite first/last two rows
    if (g == 0 || g == N - 1) {
      loadv<T, V>(R.row(i + (g == 0 ? 1 : 0)) + j, u);
      loadv<T, V>(R.row(i + (g == 0 ? 0 : -1)) + j, v);
      for (int k = 0; k < V; ++k) acc[k] = u[k] - v[k];
    } else if (g == 1 || g == N - 2) {
      loadv<T, V>(R.row(i + 1) + j, u);
      loadv<T, V>(R.row(i - 1) + j, v);
      for (int k = 0; k < V; ++k) acc[k] = (T)0.5 * (u[k] - v[k]);
    }
  } else if constexpr (OP == 7) {  // c5 rmatvec: additive
    if (g == 0) {
      loadv<T, V>(R.row(i) + j, u);
      loadv<T, V>(R.row(i + 1) + j, v);
      for (int k = 0; k < V; ++k) acc[k] -= u[k] + (T)0.5 * v[k];
    } else if (g == 1) {
      loadv<T, V>(R.row(i - 1) + j, u);
      for (int k = 0; k < V; ++k) acc[k] += u[k];
    } else if (g == 2) {
      loadv<T, V>(R.row(i - 1) + j, u);
      for (int k = 0; k < V; ++k) acc[k] += (T)0.5 * u[k];
    }
    if (g == N - 3) {
      loadv<T, V>(R.row(i + 1) + j, u);
      for (int k = 0; k < V; ++k) acc[k] -= (T)0.5 * u[k];
    } else if (g == N - 2) {
      loadv<T, V>(R.row(i + 1) + j, u);
      for (int k = 0; k < V; ++k) acc[k] -= u[k];
    } else if (g == N - 1) {
      loadv<T, V>(R.row(i - 1) + j, u);
      loadv<T, V>(R.row(i) + j, v);
      for (int k = 0; k < V; ++k) acc[k] += (T)0.5 * u[k] + v[k];
    }
  } else if constexpr (OP == 12) {  // fd2 centered matvec: overwrite
    if (g == 0) {
      loadv<T, V>(R.row(i) + j, u);
      loadv<T, V>(R.row(i + 1) + j, v);
      loadv<T, V>(R.row(i + 2) + j, w_);
      for (int k = 0; k < V; ++k) acc[k] = u[k] - (T)2 * v[k] + w_[k];
    } else if (g == N - 1) {
      loadv<T, V>(R.row(i - 2) + j, u);
      loadv<T, V>(R.row(i - 1) + j, v);
      loadv<T, V>(R.row(i) + j, w_);
      for (int k = 0; k < V; ++k) acc[k] = u[k] - (T)2 * v[k] + w_[k];
    }
  } else if constexpr (OP == 13) {  // fd2 centered rmatvec: additive
    if (g == 0) {
      loadv<T, V>(R.row(i) + j, u);
      for (int k = 0; k < V; ++k) acc[k] += u[k];
    } else if (g == 1) {
      loadv<T, V>(R.row(i - 1) + j, u);
      for (int k = 0; k < V; ++k) acc[k] -= (T)2 * u[k];
    } else if (g == 2) {
      loadv<T, V>(R.row(i - 2) + j, u);
      for (int k = 0; k < V; ++k) acc[k] += u[k];
    }
    if (g == N - 3) {
      loadv<T, V>(R.row(i + 2) + j, u);
      for (int k = 0; k < V; ++k) acc[k] += u[k];
    } else if (g == N - 2) {
      loadv<T, V>(R.row(i + 1) + j, u);
      for (int k = 0; k < V; ++k) acc[k] -= (T)2 * u[k];
    } else if (g == N - 1) {
      loadv<T, V>(R.row(i) + j, u);
      for (int k = 0; k < V; ++k) acc[k] += u[k];
    }
  }
}


#endif  // PAM_FD_DEFS_H

// CPU core for the vectorized Hungry Geese engine (host env workers).
//
// Implements the RNG-free phases of GeeseVecEnv.step() — reverse-move
// deaths, movement + food consumption, tail pop / head push, hunger
// shrink, collision deaths — as tight per-game scalar loops over the
// same struct-of-arrays numpy buffers the python engine uses (in place).
// Food REPLENISHMENT and game resets stay in python so the numpy RNG
// stream is identical in both paths: the python and native engines are
// bit-equal state machines (asserted in tests/test_vec_geese_native.py).
//
// Rules follow handyrl_amd/envs/hungry_geese.py (the single-game oracle;
// reference semantics from kaggle hungry_geese).

#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>

#include <cstdint>

namespace py = pybind11;

static constexpr int NP = 4;          // players
static constexpr int CELLS = 77;      // 7 x 11 torus
static constexpr int CAP = 77;        // body ring capacity
static constexpr int ROWS = 7, COLS = 11;
static constexpr int NFOOD = 2;
static constexpr int HUNGER_RATE = 40;

static inline int shift_cell(int cell, int a) {
    int r = cell / COLS, c = cell % COLS;
    switch (a) {
        case 0: r = (r + ROWS - 1) % ROWS; break;   // NORTH
        case 1: r = (r + 1) % ROWS; break;          // SOUTH
        case 2: c = (c + COLS - 1) % COLS; break;   // WEST
        default: c = (c + 1) % COLS; break;         // EAST
    }
    return r * COLS + c;
}

static const int OPP[4] = {1, 0, 3, 2};             // hungry_geese.OPPOSITE

void step_core(
    py::array_t<int32_t> body_a,        // (G, NP, CAP)
    py::array_t<int32_t> start_a,       // (G, NP)
    py::array_t<int32_t> length_a,      // (G, NP)
    py::array_t<bool> alive_a,          // (G, NP)
    py::array_t<int32_t> last_action_a, // (G, NP)
    py::array_t<int32_t> prev_head_a,   // (G, NP)
    py::array_t<int32_t> food_a,        // (G, NFOOD)
    py::array_t<int32_t> step_count_a,  // (G,)
    py::array_t<bool> over_a,           // (G,)
    py::array_t<uint8_t> body_grid_a,   // (G, NP, CELLS)
    py::array_t<int32_t> act_a)         // (G, NP)
{
    auto body = body_a.mutable_unchecked<3>();
    auto start = start_a.mutable_unchecked<2>();
    auto length = length_a.mutable_unchecked<2>();
    auto alive = alive_a.mutable_unchecked<2>();
    auto last_action = last_action_a.mutable_unchecked<2>();
    auto prev_head = prev_head_a.mutable_unchecked<2>();
    auto food = food_a.mutable_unchecked<2>();
    auto step_count = step_count_a.unchecked<1>();
    auto over = over_a.unchecked<1>();
    auto grid = body_grid_a.mutable_unchecked<3>();
    auto act = act_a.unchecked<2>();

    const py::ssize_t G = body_a.shape(0);
    py::gil_scoped_release release;

    for (py::ssize_t g = 0; g < G; ++g) {
        bool game_over = over(g);
        int heads[NP], new_head[NP];
        bool live[NP], ate[NP], self_crash[NP];

        // prev_head tracks ALIVE seats (finished games included, matching
        // the vectorized np.where over the full alive mask)
        for (int p = 0; p < NP; ++p) {
            int h = body(g, p, ((start(g, p) % CAP) + CAP) % CAP);
            prev_head(g, p) = alive(g, p) ? h : -1;
            heads[p] = h;
        }
        if (game_over)
            continue;

        auto kill = [&](int p) {
            alive(g, p) = false;
            length(g, p) = 0;
            for (int c = 0; c < CELLS; ++c) grid(g, p, c) = 0;
        };

        // 1) reverse-move deaths, then commit last_action for survivors
        for (int p = 0; p < NP; ++p) {
            live[p] = alive(g, p);
            int la = last_action(g, p);
            if (live[p] && la >= 0) {
                int lc = la < 0 ? 0 : (la > 3 ? 3 : la);
                if (act(g, p) == OPP[lc]) { kill(p); live[p] = false; }
            }
        }
        for (int p = 0; p < NP; ++p)
            if (live[p]) last_action(g, p) = act(g, p);

        // 2) move + food consumption (order-free: contested food implies
        // a head collision and the losers die this step anyway)
        for (int p = 0; p < NP; ++p) {
            int a = act(g, p); if (a < 0) a = 0; if (a > 3) a = 3;
            int h = heads[p]; if (h < 0) h = 0; if (h > CELLS - 1) h = CELLS - 1;
            new_head[p] = live[p] ? shift_cell(h, a) : heads[p];
            ate[p] = false;
        }
        for (int f = 0; f < NFOOD; ++f) {
            int fc = food(g, f);
            if (fc < 0) continue;
            bool any = false;
            for (int p = 0; p < NP; ++p)
                if (live[p] && new_head[p] == fc) { ate[p] = true; any = true; }
            if (any) food(g, f) = -1;
        }

        // pop tail unless fed; push new head (self-collision flagged)
        for (int p = 0; p < NP; ++p) {
            self_crash[p] = false;
            if (!live[p]) continue;
            if (!ate[p]) {
                int ti = ((start(g, p) + length(g, p) - 1) % CAP + CAP) % CAP;
                int tail = body(g, p, ti);
                length(g, p) -= 1;
                grid(g, p, tail) = 0;
            }
            int nh = new_head[p];
            self_crash[p] = grid(g, p, nh) == 1;
            int ns = ((start(g, p) - 1) % CAP + CAP) % CAP;
            start(g, p) = ns;
            body(g, p, ns) = nh;
            length(g, p) += 1;
            grid(g, p, nh) = 1;
        }

        // 3) hunger shrink every HUNGER_RATE transitions
        if ((step_count(g) + 1) % HUNGER_RATE == 0) {
            for (int p = 0; p < NP; ++p) {
                if (!live[p]) continue;
                int ti = ((start(g, p) + length(g, p) - 1) % CAP + CAP) % CAP;
                int tail = body(g, p, ti);
                length(g, p) -= 1;
                bool keep = self_crash[p] && tail == new_head[p];  // looped head
                if (!keep) grid(g, p, tail) = 0;
                if (length(g, p) <= 0) { kill(p); live[p] = false; }
            }
        }

        // 4) collision deaths: live head on a cell with >1 segment
        bool crash[NP];
        for (int p = 0; p < NP; ++p) {
            crash[p] = false;
            if (!live[p]) continue;
            int h = body(g, p, ((start(g, p) % CAP) + CAP) % CAP);
            int cnt = 0;
            for (int q = 0; q < NP; ++q) cnt += grid(g, q, h);
            crash[p] = (cnt > 1) || self_crash[p];
        }
        for (int p = 0; p < NP; ++p)
            if (crash[p]) kill(p);
    }
}

void observations_core(
    py::array_t<int32_t> body_a,        // (G, NP, CAP)
    py::array_t<int32_t> start_a,       // (G, NP)
    py::array_t<int32_t> length_a,      // (G, NP)
    py::array_t<bool> alive_a,          // (G, NP)
    py::array_t<int32_t> prev_head_a,   // (G, NP)
    py::array_t<int32_t> food_a,        // (G, NFOOD)
    py::array_t<uint8_t> body_grid_a,   // (G, NP, CELLS)
    py::array_t<uint8_t> obs_a)         // (G, 17, CELLS) canonical planes
{
    auto body = body_a.unchecked<3>();
    auto start = start_a.unchecked<2>();
    auto length = length_a.unchecked<2>();
    auto alive = alive_a.unchecked<2>();
    auto prev_head = prev_head_a.unchecked<2>();
    auto food = food_a.unchecked<2>();
    auto grid = body_grid_a.unchecked<3>();
    auto obs = obs_a.mutable_unchecked<3>();

    const py::ssize_t G = body_a.shape(0);
    py::gil_scoped_release release;

    for (py::ssize_t g = 0; g < G; ++g) {
        uint8_t* o = obs_a.mutable_data(g, 0, 0);
        // zero head/tail planes (0-7) and prev/food planes (12-16)
        for (int i = 0; i < 8 * CELLS; ++i) o[i] = 0;
        for (int i = 12 * CELLS; i < 17 * CELLS; ++i) o[i] = 0;
        for (int p = 0; p < NP; ++p) {
            if (alive(g, p)) {
                int h = body(g, p, ((start(g, p) % CAP) + CAP) % CAP);
                int ti = ((start(g, p) + length(g, p) - 1) % CAP + CAP) % CAP;
                obs(g, p, h) = 1;                       // planes 0-3: heads
                obs(g, 4 + p, body(g, p, ti)) = 1;      // planes 4-7: tails
            }
            if (prev_head(g, p) >= 0)
                obs(g, 12 + p, prev_head(g, p)) = 1;    // planes 12-15
            for (int c = 0; c < CELLS; ++c)
                obs(g, 8 + p, c) = grid(g, p, c);       // planes 8-11
        }
        for (int f = 0; f < NFOOD; ++f)
            if (food(g, f) >= 0) obs(g, 16, food(g, f)) = 1;   // plane 16
    }
}

PYBIND11_MODULE(_vec_geese_core, m) {
    m.doc() = "CPU core for the vectorized Hungry Geese engine";
    m.def("step_core", &step_core);
    m.def("observations_core", &observations_core);
}

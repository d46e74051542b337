module github.com/ashvardanian/simsimd

go 1.19
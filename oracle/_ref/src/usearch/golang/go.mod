module github.com/unum-cloud/usearch/golang

go 1.22
